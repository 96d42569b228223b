from .profile_sla import run_profile  # noqa: F401
