from byzpy_amd.engine.parameter_server.ps import ParameterServer
from byzpy_amd.engine.parameter_server.rccl import RcclParameterServer

__all__ = ["ParameterServer", "RcclParameterServer"]
