from .parser import Message, parse_line
from .watcher import Watcher
from .writer import Writer, KMSG_PATH
from .syncer import Syncer, MatchResult

__all__ = [
    "Message",
    "parse_line",
    "Watcher",
    "Writer",
    "KMSG_PATH",
    "Syncer",
    "MatchResult",
]
