CREATE TABLE le (ts TIMESTAMP TIME INDEX, name STRING PRIMARY KEY, v DOUBLE);
INSERT INTO le VALUES (1000,'alpha_1',1),(2000,'alphaX1',2),(3000,'beta%',3),(4000,'betaz',4);
SELECT name FROM le WHERE name LIKE 'alpha%' ORDER BY name;
SELECT name FROM le WHERE name LIKE 'alpha\_1' ORDER BY name;
SELECT name FROM le WHERE name LIKE 'beta\%' ORDER BY name;
SELECT name FROM le WHERE name NOT LIKE 'alpha%' ORDER BY name;
