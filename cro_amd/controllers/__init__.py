from .composableresource import ComposableResourceReconciler, ReconcileConfig  # noqa: F401
from .composabilityrequest import ComposabilityRequestReconciler  # noqa: F401
from .upstreamsyncer import UpstreamSyncer  # noqa: F401
from .setup import build_manager  # noqa: F401
