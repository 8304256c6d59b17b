from .networks import Generator, Discriminator, MappingNetwork, SynthesisNetwork  # noqa: F401
