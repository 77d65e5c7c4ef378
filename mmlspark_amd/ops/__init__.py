import torch  # noqa: F401  — loads libc10/libtorch before _hip_ops links to them
