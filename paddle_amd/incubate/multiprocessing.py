"""incubate.multiprocessing (reference: incubate/multiprocessing/ --
tensor-sharing across processes).  torch.multiprocessing provides the
same reductions over shared memory / dmabuf IPC on ROCm."""
from torch.multiprocessing import *  # noqa: F401,F403
from torch.multiprocessing import reductions  # noqa: F401
