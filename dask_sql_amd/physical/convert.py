"""RelConverter — plan-node → physical dispatch.

Mirrors the reference's physical/rel/convert.py:39-63 (dispatch keyed on
`rel.get_current_node_type()`) and utils.py Pluggable:61-91 plugin registry:
`add_plugin_class(cls, replace=True)` swaps in a converter for a node type —
the drop-in boundary of SURVEY.md §8b.
"""
from __future__ import annotations

import logging

logger = logging.getLogger(__name__)


class RelConverter:
    _plugins: dict = {}

    @classmethod
    def add_plugin_class(cls, plugin_class, replace=True):
        names = plugin_class.class_name
        if isinstance(names, str):
            names = [names]
        for name in names:
            if name in cls._plugins and not replace:
                continue
            cls._plugins[name] = plugin_class()

    @classmethod
    def convert(cls, rel, context):
        node_type = rel.get_current_node_type()
        try:
            plugin_instance = cls._plugins[node_type]
        except KeyError:
            raise NotImplementedError(
                f"No conversion for class {node_type} available (yet)"
            )
        logger.debug(
            f"Processing REL {node_type} using {plugin_instance.__class__.__name__}"
        )
        return plugin_instance.convert(rel, context=context)


class BaseRelPlugin:
    """reference physical/rel/base.py:25-124 contract."""

    class_name: str | list = None

    def convert(self, rel, context):
        raise NotImplementedError

    @staticmethod
    def assert_inputs(rel, n=1, context=None):
        # reference rel/base.py:67-86: depth-first recursion over inputs
        input_rels = rel.get_inputs()
        assert len(input_rels) == n, (len(input_rels), n)
        return [RelConverter.convert(r, context) for r in input_rels]

    @staticmethod
    def fix_column_to_row_type(cc, row_type):
        # reference rel/base.py:32-51: rename to the rel's field names
        field_names = [str(f) for f in row_type.getFieldNames()]
        cc = cc.rename(dict(zip(cc.columns, field_names)))
        return cc.limit_to(field_names)
