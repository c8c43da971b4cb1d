CREATE TABLE jm (host STRING, ts TIMESTAMP TIME INDEX, cpu DOUBLE, PRIMARY KEY (host));
CREATE TABLE jmeta (host STRING, ts TIMESTAMP TIME INDEX, team STRING, PRIMARY KEY (host));
INSERT INTO jm (host, ts, cpu) VALUES ('a', 1000, 50.0), ('b', 2000, 70.0), ('c', 3000, 90.0);
INSERT INTO jmeta (host, ts, team) VALUES ('a', 0, 'sre'), ('b', 0, 'db');
SELECT x.host, x.cpu, y.team FROM jm x JOIN jmeta y ON x.host = y.host ORDER BY x.host;
SELECT x.host, y.team FROM jm x LEFT JOIN jmeta y ON x.host = y.host ORDER BY x.host;
CREATE TABLE dv (k STRING, ts TIMESTAMP TIME INDEX, emb VECTOR(3), PRIMARY KEY (k)) WITH ('append_mode'='true');
INSERT INTO dv (k, ts, emb) VALUES ('p', 1, '[1,0,0]'), ('q', 2, '[0,1,0]'), ('r', 3, '[0.9,0.1,0]');
SELECT k, vec_l2sq_distance(emb, '[1,0,0]') AS d FROM dv ORDER BY d LIMIT 2;
