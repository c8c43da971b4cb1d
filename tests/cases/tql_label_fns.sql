CREATE TABLE tlf (ts TIMESTAMP TIME INDEX, h STRING PRIMARY KEY, v DOUBLE);
INSERT INTO tlf VALUES (30000,'web-1',5),(30000,'db-2',7);
TQL EVAL (30, 30, '30s') label_replace(tlf, 'role', '$1', 'h', '(\w+)-\d+');
TQL EVAL (30, 30, '30s') label_join(tlf, 'combo', '_', 'h', 'h');
TQL EVAL (30, 30, '30s') sum by (role) (label_replace(tlf, 'role', '$1', 'h', '(\w+)-\d+'));
