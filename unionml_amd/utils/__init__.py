from unionml_amd.utils.env import module_is_installed  # noqa: F401
