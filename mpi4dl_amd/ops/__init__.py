from .halo import TileLayout, HaloExchanger  # noqa: F401
from .spatial_conv import HaloConv2d, HaloPool2d, HaloExchangeLayer  # noqa: F401

# torchgems-compat aliases (reference src/torchgems/spatial.py)
conv_spatial = HaloConv2d
Pool = HaloPool2d
halo_exchange_layer = HaloExchangeLayer
