"""Utils: constants, names, quantities, resources, hashing, validation."""
