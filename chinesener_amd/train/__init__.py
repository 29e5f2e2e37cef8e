"""L4 training runtime: optimizers, trainer loop, metrics, checkpoints."""
