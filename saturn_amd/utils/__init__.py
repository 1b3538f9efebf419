from .ports import rendezvous_env, port_for

__all__ = ["rendezvous_env", "port_for"]
