from .pubsub import Endpoint, pub_bind, pub_connect, sub_bind, sub_connect  # noqa: F401
