from .sim import Device, Request, VLLMSim  # noqa: F401
