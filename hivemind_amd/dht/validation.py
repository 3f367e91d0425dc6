"""Pluggable DHT record validation.

Parity target: reference ``hivemind/dht/validation.py:14-122``
(``DHTRecord``, ``RecordValidatorBase`` with priority-ordered
``CompositeValidator`` and sign/strip hooks).
"""

from __future__ import annotations

import dataclasses
from typing import Iterable, List


@dataclasses.dataclass(init=True, repr=True, frozen=True)
class DHTRecord:
    key: bytes
    subkey: bytes
    value: bytes
    expiration_time: float


class DHTRecordRequestType:
    POST = "post"
    GET = "get"


class RecordValidatorBase:
    """Record validator interface: subclasses check/sign/strip stored values."""

    def validate(self, record: DHTRecord) -> bool:
        raise NotImplementedError

    def sign_value(self, record: DHTRecord) -> bytes:
        return record.value

    def strip_value(self, record: DHTRecord) -> bytes:
        return record.value

    @property
    def priority(self) -> int:
        """Validators with higher priority run earlier in a composite chain."""
        return 0

    def merge_with(self, other: "RecordValidatorBase") -> bool:
        """Try absorbing another validator of the same type; return True on success."""
        return False


class CompositeValidator(RecordValidatorBase):
    def __init__(self, validators: Iterable[RecordValidatorBase] = ()):
        self._validators: List[RecordValidatorBase] = []
        self.extend(validators)

    def extend(self, validators: Iterable[RecordValidatorBase]) -> None:
        for new_validator in validators:
            for existing in self._validators:
                if existing.merge_with(new_validator):
                    break
            else:
                self._validators.append(new_validator)
        self._validators.sort(key=lambda v: -v.priority)

    def validate(self, record: DHTRecord) -> bool:
        for i, validator in enumerate(self._validators):
            if not validator.validate(record):
                return False
            if i < len(self._validators) - 1:
                record = dataclasses.replace(record, value=validator.strip_value(record))
        return True

    def sign_value(self, record: DHTRecord) -> bytes:
        for validator in reversed(self._validators):
            record = dataclasses.replace(record, value=validator.sign_value(record))
        return record.value

    def strip_value(self, record: DHTRecord) -> bytes:
        for validator in self._validators:
            record = dataclasses.replace(record, value=validator.strip_value(record))
        return record.value

    def merge_with(self, other: RecordValidatorBase) -> bool:
        if isinstance(other, CompositeValidator):
            self.extend(other._validators)
        else:
            self.extend([other])
        return True
