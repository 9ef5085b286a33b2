from .ddp import DistributedTrainer, GradBucket  # noqa: F401
