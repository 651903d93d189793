"""modal_amd: an MI355X-native serverless function runtime.

Public API parity with the reference SDK's façade
(/root/reference/py/modal/__init__.py:12-49): App, Function/FunctionCall,
Queue, Dict, Secret, Volume, Image, Sandbox, decorators, and the dual
sync/``.aio`` calling convention — re-implemented over an in-process
scheduler that places work on the local MI355X GPUs.
"""

import sys

if sys.version_info[:2] < (3, 10):
    raise RuntimeError("modal_amd requires Python 3.10+")

__version__ = "0.1.0"

from . import billing, environments, exception, experimental, types
from ._tunnel import Tunnel, forward
from .environments import Environment
from .file_pattern_matcher import FilePatternMatcher
from .workspace import Workspace
from .app import App
from .client import Client
from .cloud_bucket_mount import CloudBucketMount
from .cls import Cls, parameter
from .config import config
from .dict import Dict
from .exception import Error
from .functions import Function, FunctionCall
from .output import enable_output
from .partial_function import (
    asgi_app,
    batched,
    concurrent,
    enter,
    exit,  # noqa: A004
    fastapi_endpoint,
    method,
    web_endpoint,
    web_server,
    wsgi_app,
)
from .image import Image
from .mount import Mount
from .network_file_system import NetworkFileSystem
from .proxy import Proxy
from .queue import Queue
from .retries import Retries
from .sandbox import ContainerProcess, FileIO, Probe, Sandbox
from .scheduler_placement import SchedulerPlacement
from .server import Server
from .snapshot import SandboxSnapshot
from .volume import FileEntry, Volume
from .runtime.execution_context import (
    current_function_call_id,
    current_input_id,
    interact,
    is_local,
)
from .schedule import Cron, Period
from .secret import Secret

__all__ = [
    "App",
    "Client",
    "CloudBucketMount",
    "Cls",
    "ContainerProcess",
    "Cron",
    "Dict",
    "Environment",
    "Error",
    "FileEntry",
    "FileIO",
    "FilePatternMatcher",
    "Function",
    "FunctionCall",
    "Image",
    "Mount",
    "NetworkFileSystem",
    "Period",
    "Probe",
    "Proxy",
    "Queue",
    "Retries",
    "Sandbox",
    "SandboxSnapshot",
    "SchedulerPlacement",
    "Secret",
    "Server",
    "billing",
    "environments",
    "Tunnel",
    "Volume",
    "Workspace",
    "experimental",
    "forward",
    "types",
    "asgi_app",
    "batched",
    "concurrent",
    "config",
    "current_function_call_id",
    "current_input_id",
    "enable_output",
    "enter",
    "exception",
    "exit",
    "fastapi_endpoint",
    "interact",
    "is_local",
    "method",
    "parameter",
    "web_endpoint",
    "web_server",
    "wsgi_app",
]
