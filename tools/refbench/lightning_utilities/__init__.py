# minimal shim of lightning_utilities for benchmarking the reference offline
from lightning_utilities.core.apply_func import apply_to_collection
from lightning_utilities.core.imports import RequirementCache, package_available
