"""Accelerator type constants for `accelerator_type=` scheduling.

Role parity: reference python/ray/util/accelerators/accelerators.py. This
build targets AMD Instinct nodes; the MI355X constant is the one that
matches real hardware here, the rest exist so configs written against the
reference import cleanly.
"""
AMD_INSTINCT_MI100 = "AMD-Instinct-MI100"
AMD_INSTINCT_MI250X = "AMD-Instinct-MI250X"
AMD_INSTINCT_MI250 = "AMD-Instinct-MI250X-MI250"
AMD_INSTINCT_MI300X = "AMD-Instinct-MI300X-OAM"
AMD_INSTINCT_MI308X = "AMD-Instinct-MI308X"
AMD_INSTINCT_MI325X = "AMD-Instinct-MI325X-OAM"
AMD_INSTINCT_MI355X = "AMD-Instinct-MI355X"
AMD_RADEON_R9_200_HD_7900 = "AMD-Radeon-R9-200-HD-7900"
AMD_RADEON_HD_7900 = "AMD-Radeon-HD-7900"
NVIDIA_TESLA_V100 = "V100"
NVIDIA_TESLA_P100 = "P100"
NVIDIA_TESLA_T4 = "T4"
NVIDIA_TESLA_P4 = "P4"
NVIDIA_TESLA_K80 = "K80"
NVIDIA_TESLA_A10G = "A10G"
NVIDIA_L4 = "L4"
NVIDIA_L40S = "L40S"
NVIDIA_A100 = "A100"
NVIDIA_H100 = "H100"
NVIDIA_H200 = "H200"
NVIDIA_B200 = "B200"
