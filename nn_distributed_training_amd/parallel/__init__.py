from .comm import Communicator, NodeLayout  # noqa: F401
