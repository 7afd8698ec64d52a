from bloombee_amd.server.server import Server  # noqa: F401
