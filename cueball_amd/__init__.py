"""cueball_amd: client-side connection pooling + DNS service discovery.

A from-scratch framework with the capabilities of node-cueball
(reference lib/index.js is the facade being mirrored here): manage a
pool of connections to a multi-node service where the nodes are listed
in DNS, with health monitoring, exponential backoff, CoDel queue-delay
shedding and rich introspection.  Built on asyncio with its own Moore
FSM runtime and DNS protocol engine.
"""

from .codel import ControlledDelay
from .connection import TcpConnection, tcp_constructor
from .connection_fsm import ClaimHandle, ConnectionSlotFSM
from .errors import (ClaimHandleMisusedError, ClaimTimeoutError,
                     ConnectionClosedError, ConnectionError_,
                     ConnectionTimeoutError, CueballError, NoBackendsError,
                     PoolFailedError, PoolStoppingError)
from .events import EventEmitter
from .fsm import FSM, FSMError
from .kang import KangServer
from .metrics import Collector, create_collector
from .pool import ConnectionPool
from .pool_monitor import monitor as pool_monitor
from .queue import Queue
from .resolver import (DNSResolver, Resolver, ResolverFSM, StaticIpResolver,
                       config_for_ip_or_domain, parse_ip_or_domain,
                       resolver_for_ip_or_domain, srv_key)
from .utils import (disable_stack_traces, enable_stack_traces,
                    stack_traces_enabled)

__version__ = "0.2.0"

__all__ = [
    "ConnectionPool",
    "ConnectionSet",
    "Resolver",
    "DNSResolver",
    "StaticIpResolver",
    "ResolverFSM",
    "resolver_for_ip_or_domain",
    "config_for_ip_or_domain",
    "parse_ip_or_domain",
    "srv_key",
    "ClaimHandle",
    "ConnectionSlotFSM",
    "ControlledDelay",
    "EventEmitter",
    "FSM",
    "FSMError",
    "Queue",
    "pool_monitor",
    "enable_stack_traces",
    "disable_stack_traces",
    "stack_traces_enabled",
    "CueballError",
    "ClaimHandleMisusedError",
    "ClaimTimeoutError",
    "NoBackendsError",
    "PoolFailedError",
    "PoolStoppingError",
    "ConnectionError_",
    "ConnectionTimeoutError",
    "ConnectionClosedError",
    "TcpConnection",
    "tcp_constructor",
    "KangServer",
    "Collector",
    "create_collector",
    "HttpAgent",
    "HttpsAgent",
]


# attach-on-demand debugging (SIGUSR2 / env toggles; the analog of the
# reference's dtrace-attach detection, lib/utils.js:59-99)
from . import debug as _debug  # noqa: E402

_debug.apply_env()
install_attach_handler = _debug.install_attach_handler
remove_attach_handler = _debug.remove_attach_handler
__all__ += ["install_attach_handler", "remove_attach_handler"]


def __getattr__(name):
    # deferred imports: the agent pulls in the HTTP client machinery
    if name in ("HttpAgent", "HttpsAgent", "PingAgent"):
        from . import agent as _agent
        return getattr(_agent, name)
    if name == "ConnectionSet":
        from .connection_set import ConnectionSet
        return ConnectionSet
    raise AttributeError("module %r has no attribute %r" % (__name__, name))
