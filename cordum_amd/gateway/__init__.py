from .app import create_app
from .auth import BasicAuthProvider, Principal, TokenBucket
