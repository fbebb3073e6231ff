from .shallow_water import ShallowWater, ModelState  # noqa: F401
