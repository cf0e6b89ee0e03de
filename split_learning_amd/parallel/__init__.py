"""Distributed split-learning runtime: transports, control plane, server, clients,
schedulers, FedAvg, clustering/selection/partition."""
