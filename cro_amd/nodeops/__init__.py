from .execs import NodeExec, LocalNodeExec, MockNodeExec, ExecError  # noqa: F401
from .kfd import GPUDevice, enumerate_gpus  # noqa: F401
from .amdgpu import AmdNodeOps, MockNodeOps, GPULoadsPresent, DriverMissing  # noqa: F401
from .cdi_spec import CDISpecWriter  # noqa: F401
