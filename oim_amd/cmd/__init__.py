"""Process entry points (reference cmd/oim-registry, cmd/oim-controller,
cmd/oim-csi-driver, cmd/oimctl)."""
