from .mamba_mixer import MambaMixer
from .selective_scan import (selective_scan, selective_scan_chunked,
                             selective_scan_ref)
