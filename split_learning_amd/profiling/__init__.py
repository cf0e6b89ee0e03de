from .layer_profiler import profile_model, network_probe, write_profiling_json  # noqa: F401
