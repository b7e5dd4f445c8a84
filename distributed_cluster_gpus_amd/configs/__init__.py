from .paper import (
    paper_scenario, single_dc_scenario, DC_GPUS_LABEL, GW_ALPHABET_LABEL,
)

__all__ = ["paper_scenario", "single_dc_scenario", "DC_GPUS_LABEL", "GW_ALPHABET_LABEL"]
