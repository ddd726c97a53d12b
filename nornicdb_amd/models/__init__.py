from .bge_m3 import BgeM3Config, BgeM3Encoder  # noqa: F401
