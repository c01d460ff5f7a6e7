from unionml_amd.utils.env import module_is_installed

if module_is_installed("bentoml"):
    from unionml_amd.services.bentoml import BentoMLService  # noqa: F401
