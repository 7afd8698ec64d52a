from bloombee_amd.models.template.generate import generate_family  # noqa: F401
