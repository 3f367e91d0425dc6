"""Schema-validated DHT records via pydantic.

Parity target: reference ``hivemind/dht/schema.py:15-185`` (``SchemaValidator``
checks stored values against a pydantic model per key prefix;
``BytesWithPublicKey`` marks subkeys that carry an ``[owner:...]`` suffix).
"""

from __future__ import annotations

import re
from typing import Any, Dict, Optional, Type

import pydantic

from ..utils.logging import get_logger
from ..utils.serializer import MSGPackSerializer
from .routing import DHTID
from .validation import DHTRecord, RecordValidatorBase

logger = get_logger(__name__)


class BytesWithPublicKey(bytes):
    """Annotation for subkeys of the form ``prefix + [owner:...]``."""

    @classmethod
    def __get_validators__(cls):
        yield cls._validate

    @classmethod
    def _validate(cls, v):
        if not isinstance(v, bytes):
            raise TypeError("bytes required")
        return v

    # pydantic v2 support
    @classmethod
    def __get_pydantic_core_schema__(cls, source_type, handler):
        from pydantic_core import core_schema

        return core_schema.no_info_plain_validator_function(cls._validate)


def conbytes(**kwargs):
    return bytes  # relaxed constraint helper for schema declarations


class SchemaValidator(RecordValidatorBase):
    """Checks that a DHT record matches a declared pydantic schema.

    Each field of the schema model corresponds to one DHT key (hashed field
    name); dictionary fields map subkey -> value type. Records for keys not in
    any schema pass through (or fail, if ``allow_extra_keys=False``).
    """

    def __init__(self, schema: Type[pydantic.BaseModel], allow_extra_keys: bool = True, prefix: Optional[str] = None):
        self._patterns_to_models: Dict[bytes, Any] = {}
        self._allow_extra_keys = allow_extra_keys
        self._add_schema(schema, prefix)

    def _add_schema(self, schema: Type[pydantic.BaseModel], prefix: Optional[str]):
        fields = schema.model_fields if hasattr(schema, "model_fields") else schema.__fields__
        for field_name, field in fields.items():
            raw_key = f"{prefix}_{field_name}" if prefix else field_name
            key_id = DHTID.generate(source=raw_key).to_bytes()
            annotation = field.annotation if hasattr(field, "annotation") else field.outer_type_
            self._patterns_to_models[key_id] = annotation

    @staticmethod
    def _is_dict_annotation(annotation) -> bool:
        return getattr(annotation, "__origin__", None) is dict

    def validate(self, record: DHTRecord) -> bool:
        model = self._patterns_to_models.get(record.key)
        if model is None:
            if self._allow_extra_keys:
                return True
            logger.debug("record key matches no schema and extra keys are disallowed")
            return False
        try:
            value = MSGPackSerializer.loads(record.value)
        except Exception:
            logger.debug("failed to deserialize record value for schema check")
            return False
        try:
            if self._is_dict_annotation(model):
                subkey_type, value_type = model.__args__
                if record.subkey:
                    try:
                        subkey = MSGPackSerializer.loads(record.subkey)
                    except Exception:
                        subkey = record.subkey
                    _check_type(subkey_type, subkey)
                _check_type(value_type, value)
            else:
                _check_type(model, value)
            return True
        except Exception as e:
            logger.debug(f"schema validation failed: {e}")
            return False

    @property
    def priority(self) -> int:
        return 5

    def merge_with(self, other: RecordValidatorBase) -> bool:
        if not isinstance(other, SchemaValidator):
            return False
        self._patterns_to_models.update(other._patterns_to_models)
        self._allow_extra_keys = self._allow_extra_keys or other._allow_extra_keys
        return True


class _CheckModelCache:
    cache: Dict[Any, Type[pydantic.BaseModel]] = {}


def _check_type(annotation, value):
    """Validate `value` against a typing annotation using a throwaway pydantic model."""
    if annotation is Any or annotation is None:
        return
    if annotation in (bytes, BytesWithPublicKey):
        if not isinstance(value, bytes):
            raise TypeError(f"expected bytes, got {type(value)}")
        return
    model = _CheckModelCache.cache.get(annotation)
    if model is None:
        model = pydantic.create_model("_Check", __config__=pydantic.ConfigDict(arbitrary_types_allowed=True), v=(annotation, ...))
        _CheckModelCache.cache[annotation] = model
    model(v=value)
