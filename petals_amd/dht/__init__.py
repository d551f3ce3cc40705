from petals_amd.dht.node import DHT, DHTNode
