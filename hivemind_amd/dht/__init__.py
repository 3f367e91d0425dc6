from .crypto import Ed25519SignatureValidator, SignatureValidator
from .dht import DHT
from .node import Blacklist, DHTNode
from .protocol import DHTProtocol
from .routing import DHTID, BinaryDHTValue, DHTKey, KBucket, RoutingTable, Subkey
from .schema import BytesWithPublicKey, SchemaValidator
from .storage import DHTLocalStorage, DictionaryDHTValue
from .traverse import simple_traverse_dht, traverse_dht
from .validation import CompositeValidator, DHTRecord, RecordValidatorBase
