"""Distributed (actor-lifecycle) callbacks (reference callback.py:14-110)."""

from typing import List, Optional, Sequence


class DistributedCallback:
    """Hooks invoked on every actor around its lifecycle events."""

    def on_init(self, actor, *args, **kwargs):
        pass

    def before_data_loading(self, actor, data, *args, **kwargs):
        pass

    def after_data_loading(self, actor, data, *args, **kwargs):
        pass

    def before_train(self, actor, *args, **kwargs):
        pass

    def after_train(self, actor, result_dict, *args, **kwargs):
        pass

    def before_predict(self, actor, *args, **kwargs):
        pass

    def after_predict(self, actor, predictions, *args, **kwargs):
        pass


class DistributedCallbackContainer:
    def __init__(self, callbacks: Optional[Sequence[DistributedCallback]]):
        self.callbacks: List[DistributedCallback] = list(callbacks or [])

    def on_init(self, actor, *args, **kwargs):
        for callback in self.callbacks:
            callback.on_init(actor, *args, **kwargs)

    def before_data_loading(self, actor, data, *args, **kwargs):
        for callback in self.callbacks:
            callback.before_data_loading(actor, data, *args, **kwargs)

    def after_data_loading(self, actor, data, *args, **kwargs):
        for callback in self.callbacks:
            callback.after_data_loading(actor, data, *args, **kwargs)

    def before_train(self, actor, *args, **kwargs):
        for callback in self.callbacks:
            callback.before_train(actor, *args, **kwargs)

    def after_train(self, actor, result_dict, *args, **kwargs):
        for callback in self.callbacks:
            callback.after_train(actor, result_dict, *args, **kwargs)

    def before_predict(self, actor, *args, **kwargs):
        for callback in self.callbacks:
            callback.before_predict(actor, *args, **kwargs)

    def after_predict(self, actor, predictions, *args, **kwargs):
        for callback in self.callbacks:
            callback.after_predict(actor, predictions, *args, **kwargs)


class EnvironmentCallback(DistributedCallback):
    """Set environment variables inside every actor process
    (reference callback.py EnvironmentCallback)."""

    def __init__(self, env_dict: dict):
        self.env_dict = dict(env_dict)

    def on_init(self, actor, *args, **kwargs):
        import os

        os.environ.update(self.env_dict)
