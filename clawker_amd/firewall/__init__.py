from .rules import EgressRulesStore, RuleKey  # noqa: F401
from .identity import IdentityAllocator  # noqa: F401
