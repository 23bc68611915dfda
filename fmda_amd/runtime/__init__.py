from .bus import MessageBus, Topic  # noqa: F401
from .streaming import FeatureRing, StreamingPredictor  # noqa: F401
from .session import MarketCalendar, MarketSession  # noqa: F401
from .producers import FeedProducers  # noqa: F401
from .assembler import StreamAssembler  # noqa: F401
