from .common import GradDivergenceProbe, Meters, build_optimizer  # noqa: F401
