from .mamba_model import MambaModel
