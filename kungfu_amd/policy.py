"""Policy framework: before/after step/epoch hooks with trained-sample
accounting.

Reference parity: srcs/python/kungfu/tensorflow/policy/{base_policy,
policy_hook}.py — policies observe training progress and drive adaptation
(batch-size scaling, cluster resize, strategy switches) from callbacks.
"""


class BasePolicy:
    def before_train(self):
        pass

    def before_epoch(self):
        pass

    def before_step(self):
        pass

    def after_step(self):
        pass

    def after_epoch(self):
        pass

    def after_train(self):
        pass


class PolicyRunner:
    """Invokes a list of policies around the training loop and accounts
    trained samples cluster-wide (reference policy_hook.py)."""

    def __init__(self, policies, batch_size):
        self.policies = list(policies)
        self.batch_size = int(batch_size)
        self.trained_samples = 0
        self.epoch = 0
        self.step = 0

    def before_train(self):
        for p in self.policies:
            p.before_train()

    def before_epoch(self):
        for p in self.policies:
            p.before_epoch()

    def before_step(self):
        for p in self.policies:
            p.before_step()

    def after_step(self):
        from kungfu_amd import size

        self.step += 1
        self.trained_samples += self.batch_size * size()
        for p in self.policies:
            p.after_step()

    def after_epoch(self):
        self.epoch += 1
        for p in self.policies:
            p.after_epoch()

    def after_train(self):
        for p in self.policies:
            p.after_train()
