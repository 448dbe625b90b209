from .seeding import set_torch_seed  # noqa: F401
