from .gps import HydraGPSConv, PerformerAttention, redraw_performer_projections
