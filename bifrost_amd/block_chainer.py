"""BlockChainer: fluent helper for chaining blocks (reference
python/bifrost/block_chainer.py surface).

    bc = bf.BlockChainer()
    bc.blocks.my_source(...)
    bc.blocks.copy('cuda')
    bc.blocks.correlate(64)
"""

import bifrost_amd.blocks as _blocks_module

__all__ = ["BlockChainer"]


class _ChainProxy(object):
    def __init__(self, chainer, module):
        self._chainer = chainer
        self._module = module

    def __getattr__(self, name):
        func = getattr(self._module, name)

        def call(*args, **kwargs):
            if self._chainer.last_block is None:
                block = func(*args, **kwargs)
            else:
                block = func(self._chainer.last_block, *args, **kwargs)
            self._chainer.last_block = block
            return block

        return call


class BlockChainer(object):
    def __init__(self):
        self.last_block = None

    @property
    def blocks(self):
        return _ChainProxy(self, _blocks_module)

    def custom(self, func):
        def call(*args, **kwargs):
            if self.last_block is None:
                block = func(*args, **kwargs)
            else:
                block = func(self.last_block, *args, **kwargs)
            self.last_block = block
            return block
        return call

    @property
    def views(self):
        import bifrost_amd.views as _views_module
        return _ChainProxy(self, _views_module)

    def print_header(self, *args, **kwargs):
        from bifrost_amd.blocks.print_header import print_header
        block = print_header(self.last_block, *args, **kwargs)
        return block
