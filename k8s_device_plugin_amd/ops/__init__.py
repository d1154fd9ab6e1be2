from .cumask import CoreMaskAllocator, cus_for_percent, hsa_cu_mask_env  # noqa: F401
