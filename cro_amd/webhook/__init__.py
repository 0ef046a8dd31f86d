from .validator import validate_composability_request, admission_validator  # noqa: F401
