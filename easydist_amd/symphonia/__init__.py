from .torch_actor import TorchDistActor, launch_actors  # noqa: F401
