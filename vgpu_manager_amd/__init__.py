"""amd-vgpu-manager: MI355X-native GPU virtualization & sharing stack
for Kubernetes (see README.md; reference parity map in SURVEY.md)."""
from .version import VERSION

__version__ = VERSION
