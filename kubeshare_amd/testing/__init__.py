"""Test fixtures the reference never had: an in-process
fake Kubernetes API server, fake GPU inventory, and CPU loopback
substrates for the native token protocol."""
